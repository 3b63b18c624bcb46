from fugue_amd.notebook.env import NotebookSetup, setup
