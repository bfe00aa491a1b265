from agentfield_amd.cli import main

main()
