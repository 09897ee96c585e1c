from .base import Action
from .actions import (CancelAction, CreateAction, DeleteAction,
                      OptimizeAction, RefreshFullAction,
                      RefreshIncrementalAction, RefreshQuickAction,
                      RestoreAction, VacuumAction, VacuumOutdatedAction)
