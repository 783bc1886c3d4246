"""Placeholder until the managed-jobs controller lands (this round)."""
def launch(task, name=None):
    raise NotImplementedError("managed jobs controller not yet wired")
def queue():
    return []
def cancel(job_ids=None, all_jobs=False):
    return 0
