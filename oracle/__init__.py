# ORACLE — TEST INFRASTRUCTURE ONLY.
# Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
# import or execute anything in this package, and only as the parity checker /
# reported CPU baseline — never as the shipped compute path. The product path
# (parseable_amd + libgpuq.so) must fail loudly when the HIP extension is
# missing; it never falls back to this code.
