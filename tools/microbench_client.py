#!/usr/bin/env python3
"""Micro-benchmark of the per-client training step on the GPU box.

Times each layer of the client hot path in isolation to locate host
overhead (bench shows ~25 ms/client for ~2 ms of GPU work):
  1. raw hipGraph replays
  2. replay + static copy-in (run_batch)
  3. dataloader iteration only
  4. full graphed epoch (run_train_epoch)
  5. full process_round
  6. eager epoch (no graphs) for comparison
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, n, sync=True):
    fn()  # warm
    if sync:
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    if sync:
        torch.cuda.synchronize()
    return (time.time() - t0) / n * 1000  # ms


def main():
    assert torch.cuda.is_available()
    from bench import build_config
    import argparse
    args = argparse.Namespace(warmup=1, steps=1, clients_per_round=10)
    config = build_config(args)
    config["model_path"] = "/tmp/mb_models"
    os.makedirs(config["model_path"], exist_ok=True)

    from msrflute_amd.core import client as client_mod
    from msrflute_amd.core.client import Client, ClientExecutor
    from msrflute_amd.models import make_model
    from msrflute_amd.models.generic_data import ArrayDataset
    from msrflute_amd.ops.arena import ParameterArena
    from tools.create_data import make_femnist_blob

    blob = make_femnist_blob(n_users=50, samples_per_user=100, seed=7)
    ds = ArrayDataset(blob, test_only=False, user_idx=-1, args={},
                      x_shape=(28, 28))
    ds.user_data = blob["user_data"]
    ds.user_data_label = blob["user_data_label"]
    client_mod.train_dataset = ds

    torch.manual_seed(1234)
    model = make_model(config["model_config"])
    server_arena = ParameterArena(model, bind_grads=True)
    ex = ClientExecutor(config, "cv_cnn_femnist", None, server_arena)

    res = {}

    # full process_round
    def full_round(i=[0]):
        c = Client([i[0] % 50], config, True)
        ex.process_round(c, 0.1, i[0])
        i[0] += 1
    res["process_round_ms"] = timeit(full_round, 30)

    # build one dataloader + graph objects directly
    from msrflute_amd.utils.dataloaders_utils import make_train_dataloader
    client_id, strcts, _, _ = Client([0], config, True).get_client_data()
    data_config = config["client_config"]["data_config"]["train"]
    dl = make_train_dataloader(data_config, None, task="cv_cnn_femnist",
                               clientx=0, data_strct=strcts[0])
    dl.to_device()

    def iterate_only():
        for b in dl:
            pass
    res["dataloader_iter_ms"] = timeit(iterate_only, 50)

    cache = ex.graph_cache
    assert cache is not None and cache.supports(), "graph path inactive!"
    batch = next(iter(dl))
    g = cache.get(batch["x"], batch["y"])

    def replays5():
        for _ in range(5):
            g.graph.replay()
    res["graph_replay5_ms"] = timeit(replays5, 50)

    def run_batch5():
        for _ in range(5):
            g.run_batch(batch["x"], batch["y"])
    res["run_batch5_ms"] = timeit(run_batch5, 50)

    def sync_loss():
        float(g.loss_acc)
    res["loss_sync_ms"] = timeit(sync_loss, 50, sync=False)

    # pure enqueue cost of a replay (no sync): is hipGraphLaunch the wall?
    res["replay_enqueue_ms"] = timeit(lambda: g.graph.replay(), 400,
                                      sync=False)
    torch.cuda.synchronize()
    from msrflute_amd.ops.graphs import epoch_graph_for
    ds_dev = dl.dataset
    eg = epoch_graph_for(cache, ds_dev.x, ds_dev.y, 20)
    if eg is not None:
        res["epoch_replay_enqueue_ms"] = timeit(
            lambda: eg.graph.replay(), 200, sync=False)
        torch.cuda.synchronize()
        order = torch.randperm(len(ds_dev.x))
        res["epoch_run_synced_ms"] = timeit(
            lambda: eg.run_epoch(ds_dev.x, ds_dev.y, order), 50)

    # graphed epoch via trainer
    from msrflute_amd.core.trainer import Trainer
    opt = ex._make_optimizer(0.1)
    tr = Trainer(model=ex.model, optimizer=opt, ss_scheduler=None,
                 train_dataloader=dl, server_replay_config=config["client_config"],
                 max_grad_norm=10.0, anneal_config=None, ignore_subtask=False,
                 arena=ex.arena)
    tr.graph_cache = cache

    def graphed_epoch():
        tr.train_desired_samples(desired_max_samples=100000)
    res["graphed_epoch_ms"] = timeit(graphed_epoch, 30)

    tr2 = Trainer(model=ex.model, optimizer=ex._make_optimizer(0.1),
                  ss_scheduler=None, train_dataloader=dl,
                  server_replay_config=config["client_config"],
                  max_grad_norm=10.0, anneal_config=None,
                  ignore_subtask=False, arena=ex.arena)
    tr2.graph_cache = None

    def eager_epoch():
        tr2.train_desired_samples(desired_max_samples=100000)
    res["eager_epoch_ms"] = timeit(eager_epoch, 30)

    # seed cost
    def reseed():
        torch.manual_seed(42)
    res["manual_seed_ms"] = timeit(reseed, 50, sync=False)

    # arena copy cost
    def copyin():
        ex.arena.copy_data_(server_arena.data)
    res["arena_copy_ms"] = timeit(copyin, 50)

    print(json.dumps(res, indent=1))


if __name__ == "__main__" and not os.environ.get("MB_THREADS"):
    main()


def thread_test():
    """Does hipGraphLaunch release the GIL? 8 epoch replays serial vs on
    4 threads."""
    import json
    from concurrent.futures import ThreadPoolExecutor
    main_res = {}
    # rebuild minimal state
    import argparse
    from bench import build_config
    from msrflute_amd.core import client as client_mod
    from msrflute_amd.core.client import Client, ClientExecutor
    from msrflute_amd.models import make_model
    from msrflute_amd.models.generic_data import ArrayDataset
    from msrflute_amd.ops.arena import ParameterArena
    from msrflute_amd.ops.graphs import epoch_graph_for
    from tools.create_data import make_femnist_blob

    args = argparse.Namespace(warmup=1, steps=1, clients_per_round=10)
    config = build_config(args)
    config["model_path"] = "/tmp/mb_models"
    os.makedirs(config["model_path"], exist_ok=True)
    blob = make_femnist_blob(n_users=8, samples_per_user=100, seed=7)
    ds = ArrayDataset(blob, test_only=False, user_idx=-1, args={},
                      x_shape=(28, 28))
    ds.user_data = blob["user_data"]
    ds.user_data_label = blob["user_data_label"]
    client_mod.train_dataset = ds

    torch.manual_seed(1)
    server_arena = ParameterArena(make_model(config["model_config"]),
                                  bind_grads=True)
    exs = [ClientExecutor(config, "cv_cnn_femnist", None, server_arena)
           for _ in range(4)]
    streams = [torch.cuda.Stream() for _ in exs]
    egs = []
    for ex in exs:
        x = torch.randn(100, 28, 28, device="cuda")
        y = torch.randint(0, 62, (100,), device="cuda")
        eg = epoch_graph_for(ex.graph_cache, x, y, 20)
        ex.graph_cache.set_lr(0.1)
        egs.append((eg, x, y))

    def one(k):
        eg, x, y = egs[k % 4]
        with torch.cuda.stream(streams[k % 4]):
            eg.run_epoch(x, y, torch.randperm(100))

    def serial8():
        for k in range(8):
            one(k)
    main_res["serial8_ms"] = timeit(serial8, 20)

    pool = ThreadPoolExecutor(max_workers=4)

    def threaded8():
        list(pool.map(one, range(8)))
    main_res["threaded8_ms"] = timeit(threaded8, 20)
    print(json.dumps(main_res))


if __name__ == "__main__" and os.environ.get("MB_THREADS"):
    thread_test()
    sys.exit(0)
