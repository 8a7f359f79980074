#!/usr/bin/env python3
"""Entry point with the reference's CLI shape: ``python main.py --feature_type ...``."""
from video_features_amd.cli import main

if __name__ == '__main__':
    main()
