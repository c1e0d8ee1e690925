"""Client identity tests (chain/identity.py).

The reference binds one ECDSA key per client
(python-sdk/bin/get_batch_accounts.sh, main.py:96); here a per-origin
HMAC key table closes the same gap: a submission claiming origin X must
carry a tag only X's key produces, so a forged origin is rejected
deterministically on every replica (VERDICT round-1 item 9).
"""
import json

import pytest

from bflc_amd.chain.client import BcosClient, CONTRACT_ADDRESS
from bflc_amd.chain.identity import (KeyTable, scores_payload,
                                     sign_with_key, update_payload)
from bflc_amd.chain.local_chain import LocalChain
from bflc_amd.chain import records
from bflc_amd.config import FLConfig


ORIGINS = [f"node_{i}" for i in range(4)]


class TestKeyTable:
    def test_sign_verify_roundtrip(self):
        kt = KeyTable(ORIGINS, seed=42)
        tag = kt.sign("update", "node_1", 7, update_payload(100, 0.25))
        assert kt.verify("update", "node_1", 7,
                         update_payload(100, 0.25), tag)

    def test_keys_are_per_origin(self):
        kt = KeyTable(ORIGINS, seed=42)
        assert kt.key("node_0") != kt.key("node_1")

    def test_wrong_key_rejected(self):
        kt = KeyTable(ORIGINS, seed=42)
        forged = sign_with_key(kt.key("node_2"), "update", "node_1", 7,
                               update_payload(100, 0.25))
        assert not kt.verify("update", "node_1", 7,
                             update_payload(100, 0.25), forged)

    def test_tampered_fields_rejected(self):
        kt = KeyTable(ORIGINS, seed=42)
        tag = kt.sign("update", "node_1", 7, update_payload(100, 0.25))
        assert not kt.verify("update", "node_1", 8,
                             update_payload(100, 0.25), tag)  # epoch
        assert not kt.verify("update", "node_1", 7,
                             update_payload(999, 0.25), tag)  # meta
        assert not kt.verify("scores", "node_1", 7,
                             update_payload(100, 0.25), tag)  # kind
        assert not kt.verify("update", "node_1", 7,
                             update_payload(100, 0.25), None)

    def test_different_seeds_different_keys(self):
        a = KeyTable(ORIGINS, seed=1)
        b = KeyTable(ORIGINS, seed=2)
        assert a.key("node_0") != b.key("node_0")

    def test_scores_payload_canonical(self):
        assert scores_payload({"b": 1.0, "a": 2.0}) == \
            scores_payload({"a": 2.0, "b": 1.0})


def make_signed_chain(tmp_path=None):
    cfg = FLConfig(client_num=4, comm_count=1, needed_update_count=2,
                   aggregate_count=1, n_features=5, n_class=2)
    keys = KeyTable([f"node_{i}" for i in range(4)], seed=cfg.seed)
    chain = LocalChain(cfg, keys=keys)
    for o in ORIGINS:
        chain.register_node(o)
    return cfg, keys, chain


def _update_json(cfg):
    dW = [[0.1] * cfg.n_class for _ in range(cfg.n_features)]
    db = [0.1] * cfg.n_class
    return records.update_record(dW, db, n_samples=10, avg_cost=0.5)


class TestSignedChain:
    def test_honest_client_accepted(self):
        cfg, keys, chain = make_signed_chain()
        c = BcosClient(chain)
        c.set_from_account_signer("node_1")  # fetches own key
        c.sendRawTransactionGetReceipt(
            CONTRACT_ADDRESS, None, "UploadLocalUpdate",
            [_update_json(cfg), 0])
        assert chain.ledger.update_count == 1

    def test_forged_origin_rejected(self):
        cfg, keys, chain = make_signed_chain()
        attacker = BcosClient(chain)
        # binds node_1's identity but holds node_3's credential
        attacker.set_from_account_signer("node_1", key=keys.key("node_3"))
        attacker.sendRawTransactionGetReceipt(
            CONTRACT_ADDRESS, None, "UploadLocalUpdate",
            [_update_json(cfg), 0])
        assert chain.ledger.update_count == 0

    def test_unsigned_rejected_when_chain_enforces(self):
        cfg, keys, chain = make_signed_chain()
        assert not chain.upload_local_update("node_1", _update_json(cfg), 0,
                                             tag=None)
        assert chain.ledger.update_count == 0

    def test_forged_scores_rejected(self):
        cfg, keys, chain = make_signed_chain()
        honest = BcosClient(chain)
        honest.set_from_account_signer("node_1")
        honest.sendRawTransactionGetReceipt(
            CONTRACT_ADDRESS, None, "UploadLocalUpdate",
            [_update_json(cfg), 0])
        scores = json.dumps({"node_1": 0.9})
        # node_0 is the committee; the attacker forges its origin
        assert not chain.upload_scores(
            "node_0", 0, scores,
            tag=sign_with_key(keys.key("node_2"), "scores", "node_0", 0,
                              scores.encode()))
        assert chain.ledger.score_count == 0

    def test_tampered_payload_rejected(self):
        cfg, keys, chain = make_signed_chain()
        up = _update_json(cfg)
        tag = sign_with_key(keys.key("node_1"), "update", "node_1", 0,
                            up.encode())
        tampered = up.replace('"n_samples": 10', '"n_samples": 9999')
        assert tampered != up
        assert not chain.upload_local_update("node_1", tampered, 0, tag=tag)
        assert chain.ledger.update_count == 0

    def test_unsigned_chain_still_works(self):
        # back-compat: a chain without a key table accepts untagged txs
        cfg = FLConfig(client_num=4, comm_count=1, needed_update_count=2,
                       aggregate_count=1)
        chain = LocalChain(cfg)
        for o in ORIGINS:
            chain.register_node(o)
        assert chain.upload_local_update("node_1", _update_json(cfg), 0)
