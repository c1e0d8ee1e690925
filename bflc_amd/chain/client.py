"""BcosClient-shaped SDK facade.

A drop-in for the surface the reference client uses
(python-sdk/main.py:13-17, 94-96, 106, 160, 198, 207, 219, 240, 245,
320): `call`, `sendRawTransactionGetReceipt`, `set_from_account_signer`,
`finish` — dispatched to a LocalChain instead of a FISCO-BCOS node over
Channel TLS. The reference main.py control flow runs unchanged in shape
against this client (see examples/run_compat_demo.py).
"""
from __future__ import annotations

from typing import Any, List, Optional, Sequence, Tuple

from bflc_amd.chain.local_chain import LocalChain

# the reference deploys the contract at this address (main.py:80)
CONTRACT_ADDRESS = "0x0000000000000000000000000000000000005006"


class BcosClient:
    """ABI-compatible local client. `contract_abi` arguments are accepted
    and ignored (dispatch is by function name, like the selector strings
    in CommitteePrecompiled.cpp:47-52)."""

    def __init__(self, chain: LocalChain) -> None:
        self._chain = chain
        self._origin: Optional[str] = None
        self._key: Optional[bytes] = None
        self._finished = False

    def _sign(self, kind: str, epoch: int, payload: str) -> Optional[bytes]:
        if self._key is None:
            return None
        from bflc_amd.chain.identity import sign_with_key
        return sign_with_key(self._key, kind, self.origin, epoch,
                             payload.encode())

    # reference main.py:96 — one chain identity per FL client. `key` is
    # the client's own HMAC credential (chain/identity.py KeyTable);
    # when the chain enforces signatures and no key is passed, the
    # client's own key is fetched from the bootstrap table (the moral
    # equivalent of reading its node_<i>.pem, get_batch_accounts.sh).
    def set_from_account_signer(self, node_id: str,
                                key: Optional[bytes] = None) -> None:
        self._origin = node_id
        if key is None and getattr(self._chain, "keys", None) is not None \
                and self._chain.keys.knows(node_id):
            key = self._chain.keys.key(node_id)
        self._key = key

    @property
    def origin(self) -> str:
        if self._origin is None:
            raise RuntimeError("no account bound: call "
                               "set_from_account_signer(node_id) first")
        return self._origin

    # ------------------------------------------------------------------
    def call(self, to_address: str, contract_abi: Any, fn_name: str,
             args: Sequence[Any] = ()) -> Tuple:
        """Read-only call (reference BcosClient.call)."""
        self._check(to_address)
        if fn_name == "QueryState":
            return self._chain.query_state(self.origin)
        if fn_name == "QueryGlobalModel":
            return self._chain.query_global_model()
        if fn_name == "QueryAllUpdates":
            return (self._chain.query_all_updates(),)
        raise ValueError(f"unknown view function {fn_name}")

    def sendRawTransactionGetReceipt(self, to_address: str,
                                     contract_abi: Any, fn_name: str,
                                     args: Sequence[Any] = ()) -> dict:
        """Signed transaction (reference BcosClient
        .sendRawTransactionGetReceipt). Returns a receipt-shaped dict."""
        self._check(to_address)
        if fn_name == "RegisterNode":
            self._chain.register_node(self.origin)
        elif fn_name == "UploadLocalUpdate":
            update, epoch = args
            self._chain.upload_local_update(
                self.origin, update, int(epoch),
                tag=self._sign("update", int(epoch), update))
        elif fn_name == "UploadScores":
            epoch, scores = args
            self._chain.upload_scores(
                self.origin, int(epoch), scores,
                tag=self._sign("scores", int(epoch), scores))
        else:
            raise ValueError(f"unknown transaction function {fn_name}")
        return {"status": "0x0", "output": "0x"}

    def finish(self) -> None:
        self._finished = True

    # ------------------------------------------------------------------
    def _check(self, to_address: str) -> None:
        if self._finished:
            raise RuntimeError("client finished")
        if to_address.lower() != CONTRACT_ADDRESS:
            raise ValueError(f"unknown contract address {to_address}")
