module github.com/agent-bom/agentbom-go

go 1.20
