"""MI355X-native model zoo.

Stands in for the reference's third-party engines (Spark MLlib classifiers,
sklearn estimators, tf.keras models — SURVEY §2.1 L4). The reference drives
engines reflectively by module path; this package is the module path the
rebuilt executor resolves, and ``translate_module_path`` maps the reference's
``tensorflow.keras...`` paths onto it so reference client scripts keep
working in shape.
"""
from typing import Optional

# tensorflow.keras module-path prefixes -> native zoo modules
_TF_TRANSLATION = {
    "tensorflow.keras.applications": "learningorchestra_amd.models.vision",
    "tensorflow.keras.models": "learningorchestra_amd.models.zoo",
    "tensorflow.keras": "learningorchestra_amd.models.zoo",
    "tensorflow": "learningorchestra_amd.models.zoo",
    "keras": "learningorchestra_amd.models.zoo",
}


def translate_module_path(module_path: str) -> Optional[str]:
    for prefix in sorted(_TF_TRANSLATION, key=len, reverse=True):
        if module_path == prefix or module_path.startswith(prefix + "."):
            return _TF_TRANSLATION[prefix]
    return None
