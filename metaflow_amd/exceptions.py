"""Exception hierarchy for metaflow_amd.

Reference parity: metaflow/exception.py (MetaflowException tree). Re-designed,
not copied: we keep only the distinctions the runtime/client actually branch on.
"""


class MFXException(Exception):
    """Base for all framework errors."""

    headline = "Flow error"

    def __init__(self, msg="", lineno=None):
        self.message = msg
        self.line_no = lineno
        super().__init__(msg)

    def __str__(self):
        prefix = "line %d: " % self.line_no if self.line_no else ""
        return "%s%s" % (prefix, self.message)


class GraphException(MFXException):
    headline = "Invalid flow graph"


class LintWarn(MFXException):
    headline = "Flow validity check failed"


class UnhandledInMergeArtifactsException(MFXException):
    headline = "Unhandled artifacts in merge"

    def __init__(self, msg, unhandled):
        super().__init__(msg)
        self.artifact_names = [k for k in unhandled]


class MissingInMergeArtifactsException(MFXException):
    headline = "Missing artifacts in merge"

    def __init__(self, msg, missing):
        super().__init__(msg)
        self.artifact_names = [k for k in missing]


class InvalidNextException(MFXException):
    headline = "Invalid self.next() transition"


class DataException(MFXException):
    headline = "Datastore error"


class DataArtifactMissingError(DataException):
    headline = "Artifact not found"


class MetadataException(MFXException):
    headline = "Metadata error"


class TaskFailedException(MFXException):
    headline = "Task failed"


class ParameterException(MFXException):
    headline = "Invalid parameter"


class NamespaceMismatchError(MFXException):
    headline = "Object not in namespace"

    def __init__(self, namespace):
        super().__init__("Object not in namespace '%s'" % namespace)


class NotFoundError(MFXException):
    headline = "Object not found"


class KernelExtensionMissing(MFXException):
    """Raised when a HIP op is requested on GPU but the native extension
    failed to load. We fail loudly instead of silently falling back to eager
    PyTorch so GPU runs always exercise the native path."""

    headline = "HIP kernel extension not loaded"
