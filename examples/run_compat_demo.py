#!/usr/bin/env python3
"""Reference-compatible BFLC demo — the python-sdk/main.py flow, TF-free.

Runs the exact reference protocol shape (reference main.py:84-358):
N client workers + 1 sponsor against the 6-function contract ABI,
each client doing  register -> poll QueryState -> local train / committee
score -> upload  with the JSON wire formats of the reference — but
against the in-process LocalChain (no FISCO-BCOS, no TLS, no solc) and
with bflc_amd's compute path instead of TensorFlow.

Differences from the reference kept deliberately:
 - workers are threads, not processes (the chain is in-process);
 - the 10-30 s poll sleeps shrink to 1-10 ms (that sleep dominated the
   reference's round wall-clock, main.py:62,231-233);
 - data is synthetic Occupancy-style tabular (no network for datasets).

Usage: python examples/run_compat_demo.py [--clients 20] [--epochs 10]
"""
import argparse
import random
import sys
import threading
import time

import numpy as np
import torch

sys.path.insert(0, __import__("os").path.dirname(
    __import__("os").path.dirname(__import__("os").path.abspath(__file__))))

from bflc_amd.chain import BcosClient, CONTRACT_ADDRESS, LocalChain, records
from bflc_amd.config import FLConfig
from bflc_amd.data import make_federated

QUERY_INTERVAL = 0.001  # reference: 10 s (main.py:62)


def local_training(client, node_index, X, Y, cfg):
    """reference main.py:103-169, numpy instead of TF1."""
    model, epoch = client.call(CONTRACT_ADDRESS, None, "QueryGlobalModel")
    model = records.deserialize(model)
    W = np.asarray(model["ser_W"], dtype=np.float32)
    b = np.asarray(model["ser_b"], dtype=np.float32)
    W0, b0 = W.copy(), b.copy()
    n = X.shape[0]
    bs = min(cfg.batch_size, n)
    lr = np.float32(cfg.learning_rate)
    avg_cost, total_batch = 0.0, max(n // bs, 1)
    for i in range(total_batch):
        xb = X[i * bs:(i + 1) * bs]
        yb = Y[i * bs:(i + 1) * bs]
        logits = xb @ W + b
        logits -= logits.max(axis=1, keepdims=True)
        e = np.exp(logits)
        p = e / e.sum(axis=1, keepdims=True)
        onehot = np.eye(cfg.n_class, dtype=np.float32)[yb]
        avg_cost += float(-np.log((p * onehot).sum(1) + 1e-12).mean()
                          ) / total_batch
        g = (p - onehot) / xb.shape[0]
        W -= lr * (xb.T @ g)
        b -= lr * g.sum(0)
    delta_W = ((W0 - W) / lr).tolist()  # main.py:153-154
    delta_b = ((b0 - b) / lr).tolist()
    update = records.update_record(delta_W, delta_b, n, avg_cost)
    client.sendRawTransactionGetReceipt(
        CONTRACT_ADDRESS, None, "UploadLocalUpdate", [update, epoch])
    return epoch


def accuracy(W, b, X, Y):
    return float(((X @ W + b).argmax(1) == Y).mean())


def local_scoring(client, X, Y, cfg):
    """reference main.py:196-229."""
    (updates,) = client.call(CONTRACT_ADDRESS, None, "QueryAllUpdates")
    if len(updates) == 0:
        return None
    updates = records.deserialize(updates)
    model, epoch = client.call(CONTRACT_ADDRESS, None, "QueryGlobalModel")
    model = records.deserialize(model)
    W0 = np.asarray(model["ser_W"], dtype=np.float32)
    b0 = np.asarray(model["ser_b"], dtype=np.float32)
    lr = np.float32(cfg.learning_rate)
    scores = {}
    for trainer_id, update in updates.items():
        up = records.parse_update(update)
        W = W0 - lr * np.asarray(up["delta_model"]["ser_W"], np.float32)
        b = b0 - lr * np.asarray(up["delta_model"]["ser_b"], np.float32)
        scores[trainer_id] = accuracy(W, b, X, Y)
    client.sendRawTransactionGetReceipt(
        CONTRACT_ADDRESS, None, "UploadScores",
        [epoch, records.serialize(scores)])
    return epoch


def run_one_node(chain, node_id, node_index, shard, cfg, max_epoch, stop):
    """reference main.py:84-277."""
    client = BcosClient(chain)
    client.set_from_account_signer(node_id)
    X = shard.x.numpy().astype(np.float32)
    Y = shard.y.numpy()
    trained_epoch = -1
    client.sendRawTransactionGetReceipt(CONTRACT_ADDRESS, None,
                                        "RegisterNode", [])
    while not stop.is_set():
        role, epoch = client.call(CONTRACT_ADDRESS, None, "QueryState")
        if epoch > max_epoch:
            break
        if epoch <= trained_epoch:
            time.sleep(random.uniform(QUERY_INTERVAL, QUERY_INTERVAL * 3))
            continue
        if role == "trainer":
            trained_epoch = local_training(client, node_index, X, Y, cfg)
        if role == "comm":
            e = local_scoring(client, X, Y, cfg)
            if e is not None:
                trained_epoch = e
        time.sleep(random.uniform(QUERY_INTERVAL, QUERY_INTERVAL * 3))
    client.finish()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--clients", type=int, default=20)
    ap.add_argument("--epochs", type=int, default=10)
    ap.add_argument("--data", default="synthetic",
                    choices=["synthetic", "occupancy"],
                    help="occupancy = the reference's real UCI CSV "
                         "(reference main.py:32-53; its published "
                         "0.9214 accuracy is on this data)")
    args = ap.parse_args()

    cfg = FLConfig() if args.clients == 20 else \
        FLConfig.for_world(args.clients)
    if args.data == "occupancy":
        from bflc_amd.data import load_occupancy
        shards, test = load_occupancy(clients=cfg.client_num)
    else:
        shards, test = make_federated(cfg)
    # per-client HMAC credentials (reference get_batch_accounts.sh):
    # the chain verifies every update/score transaction's tag
    from bflc_amd.chain.identity import KeyTable
    keys = KeyTable([f"node_{i}" for i in range(cfg.client_num)], cfg.seed)
    chain = LocalChain(cfg, log_path="bflc_chain_log.jsonl", keys=keys)

    stop = threading.Event()
    threads = []
    for i in range(cfg.client_num):
        th = threading.Thread(
            target=run_one_node,
            args=(chain, f"node_{i}", i, shards[i], cfg, args.epochs, stop),
            daemon=True)
        th.start()
        threads.append(th)

    # sponsor (reference main.py:280-340)
    sponsor = BcosClient(chain)
    sponsor.set_from_account_signer("sponsor")
    Xt = test.x.numpy().astype(np.float32)
    Yt = test.y.numpy()
    test_epoch, t0 = 0, time.time()
    while test_epoch <= args.epochs:
        model, epoch = sponsor.call(CONTRACT_ADDRESS, None,
                                    "QueryGlobalModel")
        model = records.deserialize(model)
        if epoch > test_epoch:
            W = np.asarray(model["ser_W"], np.float32)
            b = np.asarray(model["ser_b"], np.float32)
            print(f"Epoch: {test_epoch:03d}, test_acc: "
                  f"{accuracy(W, b, Xt, Yt):.4f}")  # main.py:327-328
            test_epoch = epoch
        time.sleep(QUERY_INTERVAL)
        if time.time() - t0 > 120:
            print("timeout", file=sys.stderr)
            break
    stop.set()
    for th in threads:
        th.join(timeout=5)
    print(f"done in {time.time()-t0:.2f}s "
          f"({(time.time()-t0)/max(test_epoch,1)*1e3:.0f} ms/FL-round; "
          f"the reference needs tens of seconds per round)")
    chain.close()


if __name__ == "__main__":
    main()
