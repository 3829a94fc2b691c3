#!/bin/sh
# Full Fluid113K generation driver (reference create_data.sh): 250 seeded
# scenes -> SPlisHSPlasH simulation -> 16-chunk msgpack.zst records.
# Point SPLISHSPLASH_BIN at DynamicBoundarySimulator before running
# (pass --scene-only to scene_builder.py to write scenes without it).

OUTPUT_SCENES_DIR=generate_scene
OUTPUT_DATA_DIR=generate_data
mkdir -p "$OUTPUT_SCENES_DIR" "$OUTPUT_DATA_DIR"

python scene_builder.py --output "$OUTPUT_SCENES_DIR" \
                        --seed-start 1 --num-scenes 250 \
                        --num-objects 1

python create_physics_records.py --input "$OUTPUT_SCENES_DIR" \
                                 --output "$OUTPUT_DATA_DIR"
