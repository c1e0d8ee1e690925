"""Wire-format records — key-compatible with the reference JSON schema.

The reference serializes everything as nlohmann/python JSON strings:
  Model       {"ser_W": [[...]], "ser_b": [...]}     (CommitteePrecompiled.h:24-52)
  Meta        {"n_samples": int, "avg_cost": float}  (.h:54-79)
  LocalUpdate {"delta_model": <Model json>, "meta": <Meta json>} (.h:81-107)
  scores      {trainer_id: float}                    (main.py:213-219)
Note the reference nests LocalUpdate's fields as JSON *strings* (the
`to_json_string` of each sub-struct, .h:99-104 — and main.py:155-157
builds them as objects; both appear on the wire). We accept both and
emit the object form (what python-sdk main.py:155-158 sends).
"""
from __future__ import annotations

import json
from typing import Any, Dict, List, Tuple


def serialize(data: Any) -> str:
    """reference main.py:23-26"""
    return json.dumps(data)


def deserialize(json_data: str) -> Any:
    """reference main.py:28-30"""
    return json.loads(json_data)


def model_record(ser_W: List[List[float]], ser_b: List[float]) -> str:
    return json.dumps({"ser_W": ser_W, "ser_b": ser_b})


def zero_model(n_features: int, n_class: int) -> str:
    """The on-chain initial global model (.h:30-34: zeros)."""
    return model_record([[0.0] * n_class for _ in range(n_features)],
                        [0.0] * n_class)


def parse_model(blob: str) -> Tuple[List[List[float]], List[float]]:
    d = json.loads(blob)
    return d["ser_W"], d["ser_b"]


def update_record(delta_W, delta_b, n_samples: int, avg_cost: float) -> str:
    """reference main.py:153-158"""
    return json.dumps({
        "delta_model": {"ser_W": delta_W, "ser_b": delta_b},
        "meta": {"n_samples": int(n_samples), "avg_cost": float(avg_cost)},
    })


def parse_update(blob: str) -> Dict[str, Any]:
    d = json.loads(blob)
    dm, meta = d["delta_model"], d["meta"]
    # tolerate the nested-string form of .h:99-104
    if isinstance(dm, str):
        dm = json.loads(dm)
    if isinstance(meta, str):
        meta = json.loads(meta)
    return {"delta_model": dm, "meta": meta}
