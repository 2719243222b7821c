from .docstore import DocumentStore, Collection, connect  # noqa: F401
from .metadata import Metadata  # noqa: F401
from .artifacts import ArtifactStore  # noqa: F401
from .data import Data  # noqa: F401
