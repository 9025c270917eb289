/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package commands

import (
	"github.com/spf13/cobra"

	// common imports for subcommands
	cmdgenerate "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/generate"
	cmdinit "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/init"
	cmdversion "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/version"

	// specific imports for workloads
	generateedgeplatform "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/generate/edgeplatform"
	generategateway "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/generate/gateway"
	generatemesh "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/generate/mesh"
	initedgeplatform "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/init/edgeplatform"
	initgateway "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/init/gateway"
	initmesh "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/init/mesh"
	versionedgeplatform "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/version/edgeplatform"
	versiongateway "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/version/gateway"
	versionmesh "github.com/acme/edgeplatform/cmd/edge-platform-ctl/commands/version/mesh"
	//+operator-builder:subcommands:imports
)

// EdgePlatformCtlCommand represents the base command when called without any subcommands.
type EdgePlatformCtlCommand struct {
	*cobra.Command
}

// NewEdgePlatformCtlCommand returns an instance of the EdgePlatformCtlCommand.
func NewEdgePlatformCtlCommand() *EdgePlatformCtlCommand {
	c := &EdgePlatformCtlCommand{
		Command: &cobra.Command{
			Use:   "edge-platform-ctl",
			Short: "Manage the edge platform",
			Long:  "Manage the edge platform",
		},
	}

	c.addSubCommands()

	return c
}

// Run represents the main entry point into the command
// This is called by main.main() to execute the root command.
func (c *EdgePlatformCtlCommand) Run() {
	cobra.CheckErr(c.Execute())
}

func (c *EdgePlatformCtlCommand) newInitSubCommand() {
	parentCommand := cmdinit.GetParent(cmdinit.NewBaseInitSubCommand(c.Command))
	_ = parentCommand

	// add the init subcommands
	initedgeplatform.NewEdgePlatformSubCommand(parentCommand)
	initgateway.NewEdgeGatewaySubCommand(parentCommand)
	initmesh.NewServiceMeshSubCommand(parentCommand)
	//+operator-builder:subcommands:init
}

func (c *EdgePlatformCtlCommand) newGenerateSubCommand() {
	parentCommand := cmdgenerate.GetParent(cmdgenerate.NewBaseGenerateSubCommand(c.Command))
	_ = parentCommand

	// add the generate subcommands
	generateedgeplatform.NewEdgePlatformSubCommand(parentCommand)
	generategateway.NewEdgeGatewaySubCommand(parentCommand)
	generatemesh.NewServiceMeshSubCommand(parentCommand)
	//+operator-builder:subcommands:generate
}

func (c *EdgePlatformCtlCommand) newVersionSubCommand() {
	parentCommand := cmdversion.GetParent(cmdversion.NewBaseVersionSubCommand(c.Command))
	_ = parentCommand

	// add the version subcommands
	versionedgeplatform.NewEdgePlatformSubCommand(parentCommand)
	versiongateway.NewEdgeGatewaySubCommand(parentCommand)
	versionmesh.NewServiceMeshSubCommand(parentCommand)
	//+operator-builder:subcommands:version
}

// addSubCommands adds any additional subCommands to the root command.
func (c *EdgePlatformCtlCommand) addSubCommands() {
	c.newInitSubCommand()
	c.newGenerateSubCommand()
	c.newVersionSubCommand()
}
