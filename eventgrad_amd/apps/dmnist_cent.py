"""Reference-binary-compatible entry point (dmnist-cent)."""
from ._compat import run

if __name__ == "__main__":
    raise SystemExit(run("dmnist-cent", with_trigger_args=False, with_topk=False))
