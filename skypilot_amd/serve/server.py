"""Placeholder until the serve controller lands (this round)."""
def up(task, service_name):
    raise NotImplementedError("serve controller not yet wired")
def down(service_name):
    raise NotImplementedError
def status(service_name=None):
    return []
