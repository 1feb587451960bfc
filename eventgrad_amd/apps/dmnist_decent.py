"""Reference-binary-compatible entry point (dmnist-decent)."""
from ._compat import run

if __name__ == "__main__":
    raise SystemExit(run("dmnist-decent", with_trigger_args=False, with_topk=False))
