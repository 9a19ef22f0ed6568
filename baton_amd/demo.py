"""End-to-end demo CLI — parity with /root/reference/demo.py:62-77.

Usage:
    python -m baton_amd.demo manager [--host H] [--port P]
    python -m baton_amd.demo worker  [--manager-host H] [--manager-port P] [--port P]

A manager process holds the global LinearRegressionModel; each worker
process registers, generates private synthetic data with the known true
weights (models/mlp.py), and participates in FedAvg rounds triggered via
``GET /{exp}/start_round?n_epoch=E``.
"""

from __future__ import annotations

import argparse
import logging
import random

from aiohttp import web

from baton_amd.control.manager import Manager
from baton_amd.control.worker import ExperimentWorker
from baton_amd.models.mlp import LinearRegressionModel, make_synthetic_regression
from baton_amd.utils.config import BatonConfig


class LinearTestWorker(ExperimentWorker):
    """Worker with the reference's synthetic data hook (demo.py:52-59):
    a random 5-20 batches of 32 samples per round."""

    def get_data(self):
        n = 32 * random.randint(5, 20)
        x, y = make_synthetic_regression(n, seed=random.randint(0, 2**31 - 1))
        return (x, y), n


def main() -> None:
    logging.basicConfig(level=logging.INFO)
    parser = argparse.ArgumentParser()
    parser.add_argument("role", choices=["manager", "worker"])
    parser.add_argument("--host", default="127.0.0.1")
    parser.add_argument("--port", type=int, default=8080)
    parser.add_argument("--manager-host", default="127.0.0.1")
    parser.add_argument("--manager-port", type=int, default=8080)
    args = parser.parse_args()

    config = BatonConfig()
    app = web.Application(client_max_size=1 << 30)
    model = LinearRegressionModel()

    if args.role == "manager":
        manager = Manager(app, config=config)
        manager.register_experiment(model)
    else:
        LinearTestWorker(
            app,
            model,
            manager_url=f"http://{args.manager_host}:{args.manager_port}",
            port=args.port,
            config=config,
        )
    web.run_app(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
