"""Execution / run ID formats (reference parity: utils/id_generator.go:17-29
-> exec_YYYYMMDD_HHMMSS_<8 rand>, run_...)."""
import secrets
import string
import time

_ALPHABET = string.ascii_lowercase + string.digits


def _rand(n: int = 8) -> str:
    return "".join(secrets.choice(_ALPHABET) for _ in range(n))


def _stamp() -> str:
    return time.strftime("%Y%m%d_%H%M%S")


def new_execution_id() -> str:
    return f"exec_{_stamp()}_{_rand()}"


def new_run_id() -> str:
    return f"run_{_stamp()}_{_rand()}"


def new_node_event_id() -> str:
    return f"evt_{_stamp()}_{_rand()}"
