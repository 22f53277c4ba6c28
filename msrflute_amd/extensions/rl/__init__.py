from .rl import RL, NeuralNetwork, BatchRNN

__all__ = ["RL", "NeuralNetwork", "BatchRNN"]
