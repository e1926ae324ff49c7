from .tracks import VideoStreamTrack
from .codec import select_codec, SoftwareCodec, VcnH264Codec

__all__ = ["VideoStreamTrack", "select_codec", "SoftwareCodec", "VcnH264Codec"]
