"""arroyo-amd: MI355X-native execution path for Arroyo's windowed-aggregate
operator hot path.  See DESIGN.md for the architecture and
include/arroyo_amd.h for the drop-in C ABI."""

from arroyo_amd import cabi, pipeline  # noqa: F401
