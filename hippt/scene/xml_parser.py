"""Mitsuba-like XML v1.2 scene parser (stub — implemented in milestone M5)."""
def parse_xml(path):
    raise NotImplementedError("XML scene parsing lands in a later milestone")
