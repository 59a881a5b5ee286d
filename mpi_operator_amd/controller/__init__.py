"""MPIJob v2beta1 control plane (MI355X-native rebuild of the reference Go
controller — see SURVEY.md §2.1/§3.2)."""
from .reconciler import MPIJobController, EventRecorder
from .podgroup import VolcanoCtrl, SchedulerPluginsCtrl

__all__ = ["MPIJobController", "EventRecorder", "VolcanoCtrl", "SchedulerPluginsCtrl"]
