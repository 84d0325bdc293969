#!/bin/bash
# Builds kaminpar_amd/libkaminpar_lp.so (gfx950). Run from csrc/.
set -e
cd "$(dirname "$0")"
hipcc --offload-arch=gfx950 -O3 -fPIC -std=c++17 -c lp_hip.hip -o lp_hip.o
g++ -O3 -fPIC -std=c++17 -fopenmp -Wall -c graph_gen.cpp -o graph_gen.o
g++ -O3 -fPIC -std=c++17 -fopenmp -Wall -c partition_host.cpp -o partition_host.o
hipcc -shared -fPIC lp_hip.o graph_gen.o partition_host.o -o ../libkaminpar_lp.so -lgomp -L/opt/rocm/lib -lrccl
echo "built kaminpar_amd/libkaminpar_lp.so"
