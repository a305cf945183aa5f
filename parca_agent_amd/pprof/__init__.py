from .profile import (
    DecodedProfile,
    FrameKey,
    MappingKey,
    ProfileBuilder,
    ValueType,
    decode_profile,
)

__all__ = [
    "DecodedProfile",
    "FrameKey",
    "MappingKey",
    "ProfileBuilder",
    "ValueType",
    "decode_profile",
]
