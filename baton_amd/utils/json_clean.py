"""Sanitize structures for JSON responses.

Equivalent capability to the reference's ``json_clean``
(/root/reference/utils.py:23-35): strips secrets (``key``) and tensor payloads
(``state_dict``) from dicts before they go out over HTTP, stringifies
datetimes, and recurses into containers.
"""

from __future__ import annotations

import datetime
from typing import Any

_STRIP_KEYS = frozenset({"key", "state_dict"})


def json_clean(obj: Any) -> Any:
    if isinstance(obj, dict):
        return {
            str(k): json_clean(v) for k, v in obj.items() if k not in _STRIP_KEYS
        }
    if isinstance(obj, (list, tuple, set, frozenset)):
        return [json_clean(v) for v in obj]
    if isinstance(obj, (datetime.datetime, datetime.date)):
        return obj.isoformat()
    if isinstance(obj, (str, int, float, bool)) or obj is None:
        return obj
    return str(obj)
