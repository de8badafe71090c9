from .store import Event, EventType, ObjectStore, Watch
