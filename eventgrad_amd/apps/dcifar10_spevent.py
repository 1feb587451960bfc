"""Reference-binary-compatible entry point (dcifar10-spevent)."""
from ._compat import run

if __name__ == "__main__":
    raise SystemExit(run("dcifar10-spevent", with_trigger_args=True, with_topk=True))
