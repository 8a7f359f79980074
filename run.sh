#!/usr/bin/env bash
# Example invocations (reference-parity with run.sh of the upstream repo).
set -e

# two-stream I3D with RAFT flow on 4 GPUs
python main.py --feature_type i3d --flow_type raft --device_ids 0 1 2 3 \
    --video_paths sample/v1.mp4 sample/v2.mp4 \
    --on_extraction save_numpy --output_path ./output

# CLIP features, 12 uniform frames per video, whole directory
python main.py --feature_type CLIP-ViT-B/32 --extract_method uni_12 \
    --device_ids 0 1 --video_dir ./videos --on_extraction save_pickle

# ResNet-50 per-frame features at 2 fps with big batches
python main.py --feature_type resnet50 --extraction_fps 2 --batch_size 128 \
    --device_ids 0 --video_paths sample/v1.mp4 --on_extraction save_numpy

# one long video sharded across 8 GPUs by time windows (exact)
python main.py --feature_type i3d --temporal_parallel \
    --device_ids 0 1 2 3 4 5 6 7 --video_paths long_recording.mp4 \
    --on_extraction save_numpy

# VGGish audio embeddings
python main.py --feature_type vggish --device_ids 0 \
    --video_paths sample/v1.mp4 --on_extraction print
