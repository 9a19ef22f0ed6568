from baton_amd.utils.keys import new_client_id, new_key
from baton_amd.utils.asyncio_utils import PeriodicTask, single_flight
from baton_amd.utils.progress import RunningMean
from baton_amd.utils.json_clean import json_clean

__all__ = [
    "new_client_id",
    "new_key",
    "PeriodicTask",
    "single_flight",
    "RunningMean",
    "json_clean",
]
