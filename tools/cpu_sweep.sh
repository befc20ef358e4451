#!/bin/bash
# sweep reference-CPU-baseline layouts on this host (run on the GPU box)
export MKL_THREADING_LAYER=GNU LD_LIBRARY_PATH=/opt/conda/lib
N=8192; V=512
for cfg in "4 2 2 1 64" "16 4 4 1 16" "64 8 8 1 4" "16 4 4 1 8" "8 2 2 2 32"; do
  set -- $cfg
  np=$1; px=$2; py=$3; pz=$4; omp=$5
  export OMP_NUM_THREADS=$omp
  t0=$(date +%s.%N)
  timeout 300 /opt/conda/bin/mpiexec -n $np oracle/_ref/conflux_ref $N $V $px $py $pz - /tmp/cb 1 2>/dev/null | grep _result_
  t1=$(date +%s.%N)
  echo "np=$np grid=${px}x${py}x${pz} omp=$omp wall=$(echo "$t1 $t0" | awk '{print $1-$2}')s"
done
