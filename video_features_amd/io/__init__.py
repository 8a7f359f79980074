from .sampling import (form_slices, num_samples, parse_extract_method,
                       sample_indices, timestamps_ms)
from .video import open_video
from .listing import form_list_from_user_input
from .audio import load_audio_for_video, read_wav, write_wav
from .ffmpeg import which_ffmpeg

__all__ = [
    'form_slices', 'num_samples', 'parse_extract_method', 'sample_indices',
    'timestamps_ms', 'open_video', 'form_list_from_user_input',
    'load_audio_for_video', 'read_wav', 'write_wav', 'which_ffmpeg',
]
