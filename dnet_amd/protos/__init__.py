from .wire import Frame, WireClient, WireServer, read_frame, write_frame

__all__ = ["Frame", "WireClient", "WireServer", "read_frame", "write_frame"]
