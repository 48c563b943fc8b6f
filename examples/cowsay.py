# Dependency auto-install demo: `cowsay` is not preinstalled, so the
# executor's import scan pip-installs it (from the configured wheelhouse
# / index) before running the script.
import cowsay

cowsay.cow("moo from an MI355X sandbox")
