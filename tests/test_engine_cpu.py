"""Engine CPU mode vs oracle — runs everywhere (no GPU)."""
import pytest

from layer_checks import ALL_CHECKS


@pytest.mark.parametrize("name", sorted(ALL_CHECKS))
def test_layer_cpu(name):
    ALL_CHECKS[name]("cpu")
