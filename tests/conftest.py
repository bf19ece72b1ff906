import os
import sys

import numpy as np
import pandas as pd
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X (run with -m gpu on a GPU box)")


@pytest.fixture(scope="session")
def ctx():
    from anovos_amd.shared.context import init_context

    return init_context()


@pytest.fixture(scope="session")
def income_pdf():
    """Small deterministic tabular sample mirroring the shape of the
    reference's income test dataset (numeric + categorical + nulls)."""
    rng = np.random.default_rng(42)
    n = 400
    age = rng.integers(17, 90, n).astype("float64")
    age[rng.choice(n, 20, replace=False)] = np.nan
    hours = rng.normal(40, 10, n).round(1)
    fnlwgt = rng.integers(10000, 1000000, n).astype("float64")
    wc = rng.choice(["Private", "Self-emp", "Gov", "Other"], n, p=[0.6, 0.2, 0.15, 0.05]).astype(object)
    wc[rng.choice(n, 10, replace=False)] = None
    edu = rng.choice(["HS-grad", "Bachelors", "Masters", "Doctorate", "Some-college"], n).astype(object)
    income = rng.choice(["<=50K", ">50K"], n, p=[0.75, 0.25])
    ifa = np.array([f"id_{i:05d}" for i in range(n)], dtype=object)
    return pd.DataFrame(
        {
            "ifa": ifa,
            "age": age,
            "fnlwgt": fnlwgt,
            "hours_per_week": hours,
            "workclass": wc,
            "education": edu,
            "income": income,
        }
    )


@pytest.fixture(scope="session")
def income_frame(income_pdf):
    from anovos_amd.core.frame import AnovosFrame

    return AnovosFrame.from_pandas(income_pdf)
