import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X) and the built HIP extension"
    )


@pytest.fixture(autouse=True)
def _seed_everything():
    np.random.seed(0)
    torch.manual_seed(0)


@pytest.fixture
def tiny_raw_data():
    """Three hand-built windows exercising the contract (no generator)."""
    t_register = {
        "component": "frontend", "operation": "/register", "children": [
            {"component": "user-db", "operation": "/find", "children": []},
            {"component": "user-db", "operation": "/store", "children": []},
        ],
    }
    t_compose = {
        "component": "frontend", "operation": "/compose", "children": [
            {"component": "text-svc", "operation": "/parse", "children": [
                {"component": "user-db", "operation": "/find", "children": []},
            ]},
        ],
    }
    def metrics(cpu_a, cpu_b, mem_b):
        return [
            {"component": "frontend", "resource": "cpu", "value": cpu_a},
            {"component": "user-db", "resource": "cpu", "value": cpu_b},
            {"component": "user-db", "resource": "memory", "value": mem_b},
        ]
    return [
        {"metrics": metrics(5.0, 2.0, 100.0), "traces": [t_register, t_compose]},
        {"metrics": metrics(7.5, 3.0, 110.0), "traces": [t_compose, t_compose]},
        {"metrics": metrics(1.0, 0.5, 95.0), "traces": [t_register]},
    ]
