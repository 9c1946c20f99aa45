from stoix_amd.buffers.item import ItemBuffer  # noqa: F401
from stoix_amd.buffers.trajectory import TrajectoryBuffer  # noqa: F401
from stoix_amd.buffers.per import PrioritisedBuffer  # noqa: F401
