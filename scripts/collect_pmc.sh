#!/usr/bin/env bash
# PMC counter collection recipe for round-2 kernel work (MI355X).
#
# rocprofv3 REFUSES --pmc combined with trace domains
# (-s/-r/--kernel-trace/...); collect counters in their OWN pass and
# keep kernel timing in a separate --kernel-trace --stats pass.
#
# Usage: ./scripts/collect_pmc.sh "<command>"   (run from a GPU box)
set -e
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" 2>/dev/null || cd "$(dirname "$0")/.."
mkdir -p gpurun_out/pmc
CMD=${1:-"python bench.py --steps 10 --warmup 4"}
# pass 1: timing
rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/pmc -o timing -- $CMD
# pass 2: memory + MFMA utilization counters (no trace domains!)
rocprofv3 --pmc SQ_INSTS_VALU SQ_INSTS_MFMA SQ_INSTS_LDS \
    TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum \
    --output-format csv -d gpurun_out/pmc -o pmc_mem -- $CMD
# pass 3: LDS bank conflicts
rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_BUSY_CYCLES \
    --output-format csv -d gpurun_out/pmc -o pmc_lds -- $CMD
echo "CSV outputs under gpurun_out/pmc/ — copy summaries into profiles/"
