"""``python -m video_features_amd`` — same CLI as ``python main.py``."""
from .cli import main

if __name__ == '__main__':
    main()
