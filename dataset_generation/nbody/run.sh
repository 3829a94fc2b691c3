#!/bin/sh
# Published-scale N-body generation (reference dataset_generation/nbody/run.sh):
# 5000 train trajectories of 100k isolated charged particles in 10 clusters.
# Our generator keeps the same CLI; Stick/Hinge composites via
# --n_stick/--n_hinge (rigid.py).
python -u generate_dataset.py --num-train 5000 --seed 43 \
    --n_isolated 100000 --n_workers 20 --path .. --clusters 10
