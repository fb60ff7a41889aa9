# oracle package — TEST INFRASTRUCTURE ONLY (see oracle.c header).
# Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
# import this. The product package (denormalized_amd/) must not.
