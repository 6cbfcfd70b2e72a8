#!/bin/sh
# Environment setup for an installed mlsl_amd tree (reference
# scripts/mlslvars.sh analog). Usage: source <prefix>/mlslvars.sh
MLSL_ROOT="@PREFIX@"
export MLSL_ROOT
LD_LIBRARY_PATH="${MLSL_ROOT}/lib${LD_LIBRARY_PATH:+:${LD_LIBRARY_PATH}}"
export LD_LIBRARY_PATH
PYTHONPATH="${MLSL_ROOT}/python${PYTHONPATH:+:${PYTHONPATH}}"
export PYTHONPATH
