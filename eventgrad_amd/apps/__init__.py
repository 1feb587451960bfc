"""Reference-compatible entry points (one per reference trainer binary).

Each module accepts the reference's positional CLI and maps it onto the
matching preset, e.g.

  torchrun --standalone --nproc-per-node 4 \
      -m eventgrad_amd.apps.dmnist_event 1 1 1.01

mirrors ``mpirun -np 4 ./event 1 1 1.01`` (dmnist/event/README.md:29-57).
"""
