from .settings import CHAT_MODES, FEATURES, GlobalSettings, ModelSelection, SettingsService
from .completion import (
    AutocompleteService,
    EditPredictionService,
    PredictionCache,
    build_fim_prompt,
)
from .snapshots import DiffHunk, FileSnapshot, FileSnapshotService, find_diffs
from .scm import SCMService
