"""Reference-binary-compatible entry point (dmnist-event)."""
from ._compat import run

if __name__ == "__main__":
    raise SystemExit(run("dmnist-event", with_trigger_args=True, with_topk=False))
