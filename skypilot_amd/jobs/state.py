def reconcile():
    pass
