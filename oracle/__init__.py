# ORACLE — test infrastructure only (see oracle.h).
# Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
# import this package. The product path (dgraph_amd/) must never import it.
