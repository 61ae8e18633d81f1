#!/bin/bash
# Reproduce the round's GPU evidence on an MI355X box (the commands the
# committed profiles/ artifacts came from). Run from the repo root.
set -x
python -c "import __graft_entry__ as g; g.build()"          # hipcc gfx950, in-tree .so
python -m pytest tests -m gpu -q                             # full parity suite vs oracle
python -c "import __graft_entry__ as g; g.smoke()"           # tiny reduce vs oracle
./vega_amd/host/vega_cli selftest                            # C++ host mirror goldens
python bench.py                                              # the contract line (C1, 1e9)
python bench.py --op group_count --dist zipf --steps 4 --warmup 1 --no-cpu-baseline  # C2
python bench.py --op sort  --rows 2000000000 --steps 3 --warmup 1 --no-cpu-baseline  # C3 official total size, 1 GPU
python bench.py --op join  --rows 500000000 --steps 3 --warmup 1 --no-cpu-baseline   # C4 shape
python bench.py --dtype f64 --steps 5 --warmup 2 --no-cpu-baseline                   # C1 f64 variant (deterministic)
# RCCL exchange path on hardware (1 GPU, world 1):
python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 \
  --master-port 29513 bench.py --force-dist --steps 5 --warmup 2 --no-cpu-baseline
# scatter phase breakdown (s_memtime instrumentation):
VEGA_PHASE_PROF=1 python tools/phase_prof.py --rows 200000000
# profiling (PMC in separate runs; never combined with trace domains):
export TMPDIR=/tmp
(cd /tmp && rocprofv3 --kernel-trace --stats -d $OLDPWD/gpurun_out/kt -o kt -- python $OLDPWD/bench.py --rows 200000000 --steps 2 --warmup 1 --no-cpu-baseline)
(cd /tmp && rocprofv3 --pmc FETCH_SIZE -d $OLDPWD/gpurun_out/pf -o pf -- python $OLDPWD/bench.py --rows 200000000 --steps 1 --warmup 1 --no-cpu-baseline)
(cd /tmp && rocprofv3 --pmc WRITE_SIZE -d $OLDPWD/gpurun_out/pw -o pw -- python $OLDPWD/bench.py --rows 200000000 --steps 1 --warmup 1 --no-cpu-baseline)
python tools/rocpd_summary.py gpurun_out/kt/kt_results.db
