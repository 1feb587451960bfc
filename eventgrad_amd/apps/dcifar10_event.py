"""Reference-binary-compatible entry point (dcifar10-event)."""
from ._compat import run

if __name__ == "__main__":
    raise SystemExit(run("dcifar10-event", with_trigger_args=True, with_topk=False))
