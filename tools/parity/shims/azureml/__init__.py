"""azureml namespace shim (offline) for running the reference."""
